"""`python -m pdnlp_amd` — environment / build self-check."""
import torch

import pdnlp_amd
from pdnlp_amd.ops import _try_load_ext


def main():
    print(f"pdnlp_amd {getattr(pdnlp_amd, '__version__', '0.1.0')}")
    print(f"torch {torch.__version__} | ROCm HIP {torch.version.hip}")
    ext = _try_load_ext()
    print(f"HIP extension (gfx950): {'built: ' + ext.__file__ if ext else 'NOT BUILT'}")
    if torch.cuda.is_available():
        print(f"GPUs: {torch.cuda.device_count()} x "
              f"{torch.cuda.get_device_name(0)}")
    else:
        print("GPUs: none visible (CPU/gloo test mode)")
    import torch.distributed as dist
    print(f"torch.distributed backends: nccl(RCCL)="
          f"{dist.is_nccl_available()} gloo={dist.is_gloo_available()}")


if __name__ == "__main__":
    main()
