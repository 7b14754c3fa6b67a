"""Pure-Python golden models (independent implementations used to verify the
C++ host code and the HIP kernels — SURVEY.md §5 'Unit (kernel)' tier)."""
