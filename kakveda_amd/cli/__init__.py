from kakveda_amd.cli.main import main  # noqa: F401
