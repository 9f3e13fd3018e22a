"""materialize_amd — MI355X-native incremental join/reduce engine.

A from-scratch, HIP/gfx950-native implementation of Materialize's compute
hot path (differential-dataflow join_core/reduce_core + arrangement
maintenance) behind the C ABI declared in include/mz_gpu.h, with a Python
mirror of the reference's rendering surface (render_join /
render_delta_join / render_reduce). See DESIGN.md.
"""
