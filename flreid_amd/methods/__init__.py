"""Method registry (ref:methods/__init__.py:3-14).

Each module exports Operator, Client, Server and optionally Model.
"""

import importlib

_METHOD_MODULES = {
    "baseline": "flreid_amd.methods.baseline",
    "ewc": "flreid_amd.methods.ewc",
    "mas": "flreid_amd.methods.mas",
    "icarl": "flreid_amd.methods.icarl",
    "fedavg": "flreid_amd.methods.fedavg",
    "fedprox": "flreid_amd.methods.fedprox",
    "fedcurv": "flreid_amd.methods.fedcurv",
    "fedweit": "flreid_amd.methods.fedweit",
    "fedstil": "flreid_amd.methods.fedstil",
    "fedstil-atten": "flreid_amd.methods.fedstil_atten",
}


class _LazyMethods(dict):
    """Import method modules on first access so a missing optional method
    never blocks the others."""

    def __getitem__(self, name):
        if name not in _METHOD_MODULES:
            raise KeyError(f"unknown method '{name}' "
                           f"(known: {sorted(_METHOD_MODULES)})")
        if not super().__contains__(name):
            super().__setitem__(name, importlib.import_module(_METHOD_MODULES[name]))
        return super().__getitem__(name)

    def __contains__(self, name):
        return name in _METHOD_MODULES

    def keys(self):
        return _METHOD_MODULES.keys()


methods = _LazyMethods()
