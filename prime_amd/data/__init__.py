from .loader import DataConfig, build_dataloader, SyntheticTokens, TokenFileDataset

__all__ = ["DataConfig", "build_dataloader", "SyntheticTokens", "TokenFileDataset"]
