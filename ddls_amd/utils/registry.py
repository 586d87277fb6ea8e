"""Class-path instantiation (reference ``ddls/utils.py:513-528``)."""
import importlib


def get_module_from_path(path: str):
    return importlib.import_module(path)


def get_class_from_path(path: str):
    cls_name = path.split(".")[-1]
    module_path = ".".join(path.split(".")[:-1])
    module = importlib.import_module(module_path)
    return getattr(module, cls_name)


def get_function_from_path(path: str):
    return get_class_from_path(path)
