from .local import LocalMetadataProvider


def _service_provider(flow_name, storage):
    from .service import ServiceMetadataProvider

    return ServiceMetadataProvider(flow_name, storage)


METADATA_PROVIDERS = {
    "local": LocalMetadataProvider,
    "service": _service_provider,
}

__all__ = ["LocalMetadataProvider", "METADATA_PROVIDERS"]
