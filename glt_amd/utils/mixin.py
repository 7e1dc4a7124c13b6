"""CastMixin (parity: reference python/utils/mixin.py): permissive
construction from tuples/dicts/instances."""


class CastMixin:
    @classmethod
    def cast(cls, *args, **kwargs):
        if len(args) == 1 and len(kwargs) == 0:
            x = args[0]
            if x is None:
                return None
            if isinstance(x, cls):
                return x
            if isinstance(x, (tuple, list)):
                return cls(*x)
            if isinstance(x, dict):
                return cls(**x)
        return cls(*args, **kwargs)
