"""`python -m sagecal_amd` — tool index."""
import sys

TOOLS = {
    'sagecal': 'single-node calibration (full-batch / stochastic / sim)',
    'sagecal_mpi': 'multi-band consensus calibration (torchrun ranks)',
    'buildsky': 'sky model + clusters from a FITS image (or directory)',
    'create_clusters': 'cluster file from an existing sky model',
    'restore': 'render a sky model (opt. with solutions) into FITS',
    'annotate': 'DS9 region file from sky + cluster files',
    'convert_skymodel': 'LSM <-> BBS sky-model conversion',
    'uvwriter': 'recompute MS UVW (earth or lunar frame)',
    'change_freq': 'shift an MS copy to a new centre frequency',
    'create_beam_model': 'fit element patterns into beam coefficients',
}


def main():
    if len(sys.argv) > 1 and sys.argv[1] in TOOLS:
        import importlib
        mod = importlib.import_module(f'sagecal_amd.apps.{sys.argv[1]}')
        return mod.main(sys.argv[2:])
    print("usage: python -m sagecal_amd <tool> [args]\n\ntools:")
    for k, v in TOOLS.items():
        print(f"  {k:18s} {v}")
    return 0 if len(sys.argv) == 1 else 1


if __name__ == '__main__':
    sys.exit(main())
