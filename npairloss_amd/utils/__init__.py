from .caffemodel import (
    read_caffemodel,
    write_caffemodel,
    save_caffemodel,
    load_caffemodel_into,
)

__all__ = ["read_caffemodel", "write_caffemodel", "save_caffemodel", "load_caffemodel_into"]
