"""`python -m gordo_amd` == the `gordo` CLI."""
from .cli import gordo

if __name__ == "__main__":
    gordo()
