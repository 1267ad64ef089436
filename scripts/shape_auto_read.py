"""Generate <shape> XML blocks for every OBJ in a directory.

Capability parity: reference scripts/shape_auto_read.py (XML shape-block
generation from obj dirs).

Usage: python scripts/shape_auto_read.py scene/meshes/cbox --material white
"""
import argparse
import os


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("directory")
    ap.add_argument("--material", default="white")
    ap.add_argument("--relative-to", default=None,
                    help="emit filenames relative to this dir (default: the dir itself)")
    args = ap.parse_args()
    rel = args.relative_to or os.path.dirname(args.directory.rstrip("/"))
    for fn in sorted(os.listdir(args.directory)):
        if not fn.endswith(".obj"):
            continue
        path = os.path.relpath(os.path.join(args.directory, fn), rel)
        print(f'    <shape type="obj">\n'
              f'        <string name="filename" value="{path}"/>\n'
              f'        <ref type="material" id="{args.material}"/>\n'
              f'    </shape>')


if __name__ == "__main__":
    main()
