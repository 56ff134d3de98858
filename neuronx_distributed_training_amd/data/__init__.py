from .datamodule import (  # noqa: F401
    BaseDataModule,
    HFDataModule,
    SyntheticDataModule,
    build_datamodule,
)
