"""CDNA4 HIP kernel layer (MI355X/gfx950).

`kernels` — GPU launch wrappers (fail loudly without the extension);
`reference` — plain-PyTorch fp32 oracle used by parity tests;
`_build` — hipcc build driver producing the in-tree libinfomesh_hip.so.
"""
from . import _ext

def extension_available() -> bool:
    return _ext.available()
