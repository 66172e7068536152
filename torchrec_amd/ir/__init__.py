"""IR export (reference: torchrec/ir/__init__.py)."""

from torchrec_amd.ir.serializer import (  # noqa: F401
    decapsulate_ir_modules,
    encapsulate_ir_modules,
)
