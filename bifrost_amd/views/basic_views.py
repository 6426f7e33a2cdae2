"""Basic header-transform views (reference views/basic_views.py
semantics): each returns a shallow block copy whose output rings apply
a header transform on the fly (pipeline.block_view)."""

import numpy as np

from bifrost_amd.DataType import DataType
from bifrost_amd.pipeline import block_view
from bifrost_amd.units import convert_units

__all__ = ["custom", "rename_axis", "reinterpret_axis", "reverse_scale",
           "add_axis", "delete_axis", "astype", "split_axis",
           "merge_axes"]


def custom(block, hdr_transform):
    """Alias for `bifrost_amd.pipeline.block_view`."""
    return block_view(block, hdr_transform)


def rename_axis(block, old, new):
    def header_transform(hdr, old=old, new=new):
        axis = hdr["_tensor"]["labels"].index(old)
        hdr["_tensor"]["labels"][axis] = new
        return hdr
    return block_view(block, header_transform)


def reinterpret_axis(block, axis, label, scale=None, units=None):
    """Manually reinterpret the label/scale/units of an axis."""
    def header_transform(hdr, axis=axis, label=label, scale=scale,
                         units=units):
        tensor = hdr["_tensor"]
        if isinstance(axis, str):
            axis = tensor["labels"].index(axis)
        if label is not None:
            tensor["labels"][axis] = label
        if scale is not None:
            # copy: the header may be transformed repeatedly and later
            # transforms mutate scale lists in place
            tensor["scales"][axis] = list(scale)
        if units is not None:
            tensor["units"][axis] = units
        return hdr
    return block_view(block, header_transform)


def reverse_scale(block, axis):
    """Manually reverse the scale step of an axis."""
    def header_transform(hdr, axis=axis):
        tensor = hdr["_tensor"]
        if isinstance(axis, str):
            axis = tensor["labels"].index(axis)
            tensor["scales"][axis][1] *= -1
        return hdr
    return block_view(block, header_transform)


def add_axis(block, axis, label=None, scale=None, units=None):
    """Insert a length-1 dimension at `axis` (after the named axis when
    a string is given)."""
    def header_transform(hdr, axis=axis, label=label, scale=scale,
                         units=units):
        tensor = hdr["_tensor"]
        if isinstance(axis, str):
            axis = tensor["labels"].index(axis) + 1
        if axis < 0:
            axis += len(tensor["shape"]) + 1
        tensor["shape"].insert(axis, 1)
        if "labels" in tensor:
            tensor["labels"].insert(axis, label)
        if "scales" in tensor:
            tensor["scales"].insert(axis,
                                    list(scale) if scale is not None
                                    else None)
        if "units" in tensor:
            tensor["units"].insert(axis, units)
        return hdr
    return block_view(block, header_transform)


def delete_axis(block, axis):
    """Remove a length-1 dimension at `axis`."""
    def header_transform(hdr, axis=axis):
        tensor = hdr["_tensor"]
        specified = axis
        if isinstance(axis, str):
            axis = tensor["labels"].index(axis)
        if axis < 0:
            axis += len(tensor["shape"]) + 1
        if tensor["shape"][axis] != 1:
            raise ValueError("Cannot delete non-unitary axis %r with "
                             "shape %s" % (specified,
                                           tensor["shape"][axis]))
        del tensor["shape"][axis]
        for key in ("labels", "scales", "units"):
            if key in tensor:
                del tensor[key][axis]
        return hdr
    return block_view(block, header_transform)


def astype(block, dtype):
    """Reinterpret the data type (adjusting the fastest dim size)."""
    def header_transform(hdr, new_dtype=dtype):
        tensor = hdr["_tensor"]
        old_itemsize = DataType(tensor["dtype"]).itemsize
        new_itemsize = DataType(new_dtype).itemsize
        old_axissize = old_itemsize * tensor["shape"][-1]
        if old_axissize % new_itemsize:
            raise ValueError("New type not compatible with data shape")
        tensor["shape"][-1] = old_axissize // new_itemsize
        tensor["dtype"] = str(DataType(new_dtype))
        return hdr
    return block_view(block, header_transform)


def split_axis(block, axis, n, label=None):
    """Split `axis` into (axis, n) pairs of dims."""
    def header_transform(hdr, axis=axis, n=n, label=label):
        tensor = hdr["_tensor"]
        if isinstance(axis, str):
            axis = tensor["labels"].index(axis)
        shape = tensor["shape"]
        if shape[axis] == -1:  # frame axis
            hdr["gulp_nframe"] = (hdr["gulp_nframe"] - 1) // n + 1
        else:
            if shape[axis] % n:
                raise ValueError("Split does not evenly divide axis "
                                 "(%s // %s)" % (shape[axis], n))
            shape[axis] //= n
        shape.insert(axis + 1, n)
        if "units" in tensor:
            tensor["units"].insert(axis + 1, tensor["units"][axis])
        if "labels" in tensor:
            if label is None:
                label = tensor["labels"][axis] + "_split"
            tensor["labels"].insert(axis + 1, label)
        if "scales" in tensor:
            tensor["scales"].insert(axis + 1,
                                    [0, tensor["scales"][axis][1]])
            tensor["scales"][axis][1] *= n
        return hdr
    return block_view(block, header_transform)


def merge_axes(block, axis1, axis2, label=None):
    """Merge two adjacent dims into one."""
    def header_transform(hdr, axis1=axis1, axis2=axis2, label=label):
        tensor = hdr["_tensor"]
        if isinstance(axis1, str):
            axis1 = tensor["labels"].index(axis1)
        if isinstance(axis2, str):
            axis2 = tensor["labels"].index(axis2)
        axis1, axis2 = sorted([axis1, axis2])
        if axis2 != axis1 + 1:
            raise ValueError("Merge axes must be adjacent")
        n = tensor["shape"][axis2]
        if n == -1:
            raise ValueError("Second merge axis cannot be frame axis")
        elif tensor["shape"][axis1] == -1:
            hdr["gulp_nframe"] *= n
        else:
            tensor["shape"][axis1] *= n
        del tensor["shape"][axis2]
        if "scales" in tensor and "units" in tensor:
            scale1 = tensor["scales"][axis1][1]
            scale2 = tensor["scales"][axis2][1]
            scale2 = convert_units(scale2, tensor["units"][axis2],
                                   tensor["units"][axis1])
            if not np.isclose(scale1, n * scale2):
                raise ValueError("Scales of merge axes do not line up: "
                                 "%s != %s" % (scale1, n * scale2))
            tensor["scales"][axis1][1] = scale2
            del tensor["scales"][axis2]
            del tensor["units"][axis2]
        if "labels" in tensor:
            if label is not None:
                tensor["labels"][axis1] = label
            del tensor["labels"][axis2]
        return hdr
    return block_view(block, header_transform)
