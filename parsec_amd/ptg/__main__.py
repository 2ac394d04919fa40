"""`python -m parsec_amd.ptg <file.jdf>` — the parsec_ptgpp driver.

Counterpart of the reference's ptg-compiler/main.c CLI: compiles a .jdf
into a loadable gfx950 module (cached by content hash) and prints the
paths. Options:
  --emit-cpp   print the generated C++ to stdout instead of compiling
  -v           verbose
"""
import argparse
import sys


def main():
    ap = argparse.ArgumentParser(prog="parsec_ptgpp")
    ap.add_argument("jdf", help=".jdf source file")
    ap.add_argument("--emit-cpp", action="store_true",
                    help="print generated C++ and exit")
    ap.add_argument("-v", "--verbose", action="store_true")
    args = ap.parse_args()

    from parsec_amd.ptg import compile_jdf, generate_cpp, parse_jdf
    if args.emit_cpp:
        import os
        import re
        with open(args.jdf) as f:
            text = f.read()
        name = re.sub(r"\W", "_",
                      os.path.splitext(os.path.basename(args.jdf))[0])
        sys.stdout.write(generate_cpp(parse_jdf(text), name))
        return
    mod = compile_jdf(args.jdf, verbose=args.verbose)
    print(mod.so_path)


if __name__ == "__main__":
    main()
