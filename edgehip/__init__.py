"""edgehip — MI355X-native FUSE object-storage data path.

The product is native (C++/HIP): the `edgehip` CLI mounts the filesystem and
libedgehip.so carries the data path. This Python package is the test/bench
harness only (ctypes bindings + golden models + mount control); nothing in the
data path runs in Python.
"""

from .lib import EhLib, get_lib  # noqa: F401
from .mount import Mount  # noqa: F401
