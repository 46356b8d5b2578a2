from ._operation import all_to_all_uneven

__all__ = ["all_to_all_uneven"]
