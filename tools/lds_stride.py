import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch
from agentfield_amd.ops import _lib

src = (torch.arange(256, dtype=torch.uint8, device="cuda") + 1)
for size in (1, 2, 4):
    out = torch.zeros(1024, dtype=torch.uint8, device="cuda")
    _lib.check(_lib.lib().af_lds_stride_probe(
        _lib.ptr(out), _lib.ptr(src), size, _lib.cur_stream()), "probe")
    torch.cuda.synchronize()
    o = out.cpu()
    nz = (o != 0xEE).nonzero().flatten()
    print(f"size={size}: wrote {len(nz)} bytes; first 16 idx/val:",
          [(int(i), int(o[i])) for i in nz[:16]])
