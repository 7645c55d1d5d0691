"""``python -m clearml_serving_amd`` == the clearml-serving-amd CLI."""

from .cli import main

if __name__ == "__main__":
    main()
