from kungfu_amd.launcher.run import main as kungfu_run_main  # noqa
