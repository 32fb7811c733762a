from .mnist_data import (DataSet, Datasets, SyntheticDataSet, load_mnist,  # noqa: F401
                         read_data_sets)
