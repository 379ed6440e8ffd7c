from .training import RegressionDataset, RegressionModel
