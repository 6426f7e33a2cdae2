"""Header-transform views (reference python/bifrost/views/basic_views.py
surface): wrap a block so downstream blocks see a transformed sequence
header — no data movement."""

from bifrost_amd.views.basic_views import (astype, split_axis,  # noqa: F401
                                           merge_axes, rename_axis,
                                           delete_axis, add_axis, custom,
                                           reinterpret_axis, reverse_scale)
