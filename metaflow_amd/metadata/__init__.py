from .local import LocalMetadataProvider, METADATA_PROVIDERS

__all__ = ["LocalMetadataProvider", "METADATA_PROVIDERS"]
