"""`python -m kungfu_amd.run` == kungfu-run."""
from kungfu_amd.launcher.run import main

if __name__ == "__main__":
    main()
