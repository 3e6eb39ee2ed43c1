"""Batch-element containers (reference: libai/data/structures.py:32-196).

``DistTensorData`` keeps the reference's metadata contract — which pipeline
stage a field belongs to (placement_idx: 0 = first, -1 = last) — but in the
explicit-mesh world it is just a tagged tensor: every rank loads the batch
and the pipeline engine reads only the fields its stage needs.
"""

import torch

__all__ = ["DistTensorData", "Instance"]


class DistTensorData:
    def __init__(self, tensor, sbp_list=None, placement_idx=0):
        self.tensor = tensor
        self.sbp_list = sbp_list or ["split_batch"]
        self.placement_idx = placement_idx

    def to(self, *args, **kwargs):
        self.tensor = self.tensor.to(*args, **kwargs)
        return self

    @property
    def shape(self):
        return self.tensor.shape

    def __repr__(self):
        return f"DistTensorData(shape={tuple(self.tensor.shape)}, placement_idx={self.placement_idx})"

    @staticmethod
    def stack(items):
        assert all(isinstance(x, DistTensorData) for x in items)
        t = torch.stack([x.tensor for x in items])
        return DistTensorData(t, items[0].sbp_list, items[0].placement_idx)


class Instance:
    """Named-field sample container with batch ``stack`` (reference:
    structures.py:108-196)."""

    def __init__(self, **kwargs):
        self._fields = {}
        for k, v in kwargs.items():
            self.set(k, v)

    def set(self, name, value):
        self._fields[name] = value

    def get(self, name):
        return self._fields[name]

    def has(self, name):
        return name in self._fields

    def remove(self, name):
        del self._fields[name]

    def get_fields(self):
        return self._fields

    def __setattr__(self, name, val):
        if name.startswith("_"):
            super().__setattr__(name, val)
        else:
            self.set(name, val)

    def __getattr__(self, name):
        if name.startswith("_"):
            raise AttributeError(name)
        try:
            return self._fields[name]
        except KeyError:
            raise AttributeError(f"Instance has no field {name!r}") from None

    def __contains__(self, name):
        return name in self._fields

    def __len__(self):
        return len(self._fields)

    def __repr__(self):
        return f"Instance(fields={list(self._fields.keys())})"

    @staticmethod
    def stack(instances):
        """Stack a list of Instances into one batched Instance."""
        assert len(instances) > 0
        batched = Instance()
        for key in instances[0].get_fields():
            values = [inst.get(key) for inst in instances]
            v0 = values[0]
            if isinstance(v0, DistTensorData):
                batched.set(key, DistTensorData.stack(values))
            elif torch.is_tensor(v0):
                batched.set(key, torch.stack(values))
            else:
                batched.set(key, values)
        return batched

    def to_dict(self):
        """Tensors-by-name view for model(**batch) calls."""
        out = {}
        for k, v in self._fields.items():
            out[k] = v.tensor if isinstance(v, DistTensorData) else v
        return out
