"""`python -m hippt scene.xml` == the offline render CLI (reference `pt`)."""
from .cli import main

if __name__ == "__main__":
    main()
