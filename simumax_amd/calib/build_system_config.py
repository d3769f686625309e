"""One-click MI355X system-config builder.

Parity target: tools/b200/build_current_machine_system_config.py: run the
compute sweeps (GEMM / grouped / SDP / bandwidth) on this GPU, fold into
configs/system/mi355x.json, and — when launched under torchrun with >1
rank — the RCCL collective sweeps too.

Usage (single GPU):   python -m simumax_amd.calib.build_system_config
Usage (8-GPU node):   python -m torch.distributed.run --standalone \
                        --nproc-per-node 8 -m simumax_amd.calib.build_system_config
"""

import os
import sys


def main():
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if world > 1:
        from . import rccl_sweep

        rccl_sweep.main()
        if rank != 0:
            return
    if rank == 0:
        from . import sweeps

        sys.argv = ["sweeps", "all"]
        sweeps.main()
        from . import merge

        merge.main()
        if world > 1:
            from . import merge_rccl

            merge_rccl.main()


if __name__ == "__main__":
    main()
