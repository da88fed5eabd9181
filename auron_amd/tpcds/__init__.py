from .schema import SCHEMAS, table_schema  # noqa: F401
from .datagen import generate_table, write_dataset, dataset_paths  # noqa: F401
from .queries import QUERIES  # noqa: F401
