from .ledger import InvocationLedger, InvocationRecord  # noqa: F401
