from rocps.checkpoint.bundle import (  # noqa: F401
    write_bundle, read_bundle, list_bundle, BundleWriter, BundleReader,
    TF_DTYPE, TORCH_DTYPE)
