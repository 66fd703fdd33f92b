"""Diagnostics: ``python -m torchstore_amd`` prints the environment the
transport auto-selection will see on this machine."""

import os
import socket

import torch


def main() -> None:
    import torchstore_amd
    from torchstore_amd.ops import gpu
    from torchstore_amd.transport.base import _env_on

    print(f"torchstore_amd {torchstore_amd.__version__}")
    print(f"  host: {os.environ.get('HOSTNAME') or socket.gethostname()}")
    print(f"  torch {torch.__version__} (hip {torch.version.hip})")
    print(f"  gpus visible: {torch.cuda.device_count() if torch.cuda.is_available() else 0}")
    try:
        ext = gpu.ext()
        print(f"  _hipstore extension: loaded ({ext.__file__})")
        print(f"  hip devices (ext): {ext.device_count()}")
    except Exception as exc:  # noqa: BLE001
        print(f"  _hipstore extension: NOT available ({exc})")
    gates = {
        "HIP_IPC": _env_on("TORCHSTORE_AMD_IPC_ENABLED"),
        "SHARED_MEMORY": _env_on("TORCHSTORE_AMD_SHM_ENABLED"),
        "RCCL": _env_on("TORCHSTORE_AMD_RCCL_ENABLED"),
        "GLOO": _env_on("TORCHSTORE_AMD_GLOO_ENABLED"),
    }
    print("  transport gates:", ", ".join(
        f"{k}={'on' if v else 'OFF'}" for k, v in gates.items()
    ))
    chunk_mb = os.environ.get("TORCHSTORE_AMD_IPC_CHUNK_MB", "512")
    print(f"  ipc staging chunk: {chunk_mb} MB; "
          f"sd pipeline: {os.environ.get('TORCHSTORE_AMD_SD_PIPELINE', '8')}; "
          f"mutable shm: {os.environ.get('TORCHSTORE_AMD_MUTABLE_SHM', '0')}")
    if torch.cuda.is_available():
        free, cap = torch.cuda.mem_get_info()
        print(f"  HBM: {free / 1e9:.0f} GB free / {cap / 1e9:.0f} GB")


if __name__ == "__main__":
    main()
