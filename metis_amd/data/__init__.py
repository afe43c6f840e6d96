from metis_amd.data.dataset import TokenDataset, TokenLoader  # noqa: F401
