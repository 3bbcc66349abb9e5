"""Restricted (de)serialization for control-plane RPC.

The reference pickles message dataclasses and guards unpickling with an
allow-list (ref: dlrover/python/util/dlrover_pickle.py,
dlrover/python/common/comm.py:105). We do the same: only classes from
modules on the allow-list may be instantiated from the wire.
"""

import io
import pickle

_ALLOWED_MODULE_PREFIXES = (
    "dlrover_amd.common.comm",
    "dlrover_amd.common.node",
    "dlrover_amd.diagnosis",
    "collections",
    "datetime",
)

# builtins must be NAME-allowlisted, not module-allowlisted: a blanket
# "builtins" entry would let a REDUCE opcode resolve builtins.eval/exec/
# getattr — remote code execution from the control-plane wire
_SAFE_BUILTINS = frozenset(
    {
        "set",
        "frozenset",
        "complex",
        "bytearray",
        "bytes",
        "list",
        "tuple",
        "dict",
        "int",
        "float",
        "bool",
        "str",
        "slice",
        "range",
    }
)


class _RestrictedUnpickler(pickle.Unpickler):
    def find_class(self, module, name):
        if module == "builtins":
            if name in _SAFE_BUILTINS:
                return super().find_class(module, name)
            raise pickle.UnpicklingError(
                f"dlrover_amd RPC refuses builtins.{name}"
            )
        if any(module == p or module.startswith(p + ".") for p in _ALLOWED_MODULE_PREFIXES):
            return super().find_class(module, name)
        raise pickle.UnpicklingError(
            f"dlrover_amd RPC refuses to unpickle {module}.{name}: "
            "module not on the control-plane allow-list"
        )


def dumps(obj) -> bytes:
    return pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)


def loads(data: bytes):
    return _RestrictedUnpickler(io.BytesIO(data)).load()
