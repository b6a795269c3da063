"""FunctionTransformer helpers (spec: gordo/machine/model/transformer_funcs/general.py).

Example config use:
    sklearn.preprocessing.FunctionTransformer:
        func: gordo_amd.machine.model.transformer_funcs.general.multiply_by
        kw_args:
            factor: 2
"""


def multiply_by(X, factor):
    """
    >>> multiply_by(2, 3)
    6
    """
    return X * factor
