"""Empirically map ds_read_tr16_b64: which LDS element reaches which
lane/slot, for several address patterns."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from torch.utils import cpp_extension

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
os.makedirs("/tmp/tr_probe_build", exist_ok=True)
mod = cpp_extension.load(
    name="tr_probe_ext",
    sources=[os.path.join(os.path.dirname(__file__), "..", "ray_amd",
                          "csrc", "hip", "tr_probe.hip")],
    build_directory="/tmp/tr_probe_build",
    verbose=False,
)

def run(label, addr_fn):
    addrs = torch.tensor([addr_fn(l) for l in range(64)],
                         dtype=torch.int32, device="cuda")
    out = mod.tr_probe(addrs).cpu().view(64, 4)
    print(f"== {label}")
    for l in range(0, 64, 1):
        print(f"lane {l:2d} addr {int(addrs[l]):4d} -> "
              + " ".join(f"{int(x):4d}" for x in out[l]))

run("m156: (l&15) + (l>>4)*64", lambda l: (l & 15) + (l >> 4) * 64)
run("linear: 4*l", lambda l: 4 * l)
