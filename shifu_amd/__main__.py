"""`python -m shifu_amd` — alias for the training CLI (shifu_amd.run)."""
from shifu_amd.run import main

if __name__ == "__main__":
    raise SystemExit(main())
