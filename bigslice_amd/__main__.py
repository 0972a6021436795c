"""`python -m bigslice_amd`: environment and build info."""


import torch

import bigslice_amd as bs
from bigslice_amd import kernels


def main():
    print(f"bigslice_amd {bs.__version__}")
    print(f"torch {torch.__version__} (hip {torch.version.hip})")
    print(f"HIP extension loaded: {kernels.have_extension()}")
    print(f"GPU available: {torch.cuda.is_available()}")
    if torch.cuda.is_available():
        print(f"device: {torch.cuda.get_device_name(0)}")
    print(f"registered funcs: {len(bs.runtime.session._funcs)}")


if __name__ == "__main__":
    main()
