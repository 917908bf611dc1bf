from .pyfunc import PythonModel, PyFuncModel, log_model, load_model, predict_udf, predict_table

__all__ = ["PythonModel", "PyFuncModel", "log_model", "load_model", "predict_udf", "predict_table"]
