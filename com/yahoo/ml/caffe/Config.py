"""Config facade (reference python Config.py: attribute get/set proxied
to the Scala Config via `name_$eq`; here it wraps the native Config
directly with the same constructor shape `Config(sc, args)`)."""

from caffeonspark_amd.api.config import Config as _NativeConfig


class Config:
    def __init__(self, sc=None, args=None):
        self.sc = sc
        self._native = _NativeConfig(list(args or []))

    def __getattr__(self, name):
        if name.startswith("_") or name == "sc":
            raise AttributeError(name)
        return getattr(self._native, name)

    def __setattr__(self, name, value):
        if name in ("sc", "_native"):
            object.__setattr__(self, name, value)
        else:
            setattr(self._native, name, value)

    @property
    def native(self):
        return self._native
