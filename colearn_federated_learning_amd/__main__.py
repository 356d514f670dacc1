"""``python -m colearn_federated_learning_amd`` == the cli entry point."""

from .cli import main

main()
