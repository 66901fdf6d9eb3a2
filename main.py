"""Drop-in root entrypoint matching the reference layout
(/root/reference/main.py): `python main.py [flags]` from the repo root
trains exactly as `python -m factorvae_amd.main [flags]` does.
"""
from factorvae_amd.main import build_argparser, main

if __name__ == "__main__":
    main()
