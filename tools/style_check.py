"""Style gate (reference tools/style_check.py:22-26 ran pycodestyle over
the core, excluding examples).  pycodestyle is not available offline, so
this enforces the load-bearing subset directly: line length <= 79 for
core files (<= 100 for tests/scripts), no trailing whitespace, no tabs,
and every core module compiles."""

import os
import py_compile
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CORE = os.path.join(REPO, 'parallax_amd')


def check_file(path, max_len):
    errors = []
    with open(path) as f:
        for i, line in enumerate(f, 1):
            raw = line.rstrip('\n')
            if len(raw) > max_len:
                errors.append('%s:%d: line too long (%d > %d)'
                              % (path, i, len(raw), max_len))
            if raw != raw.rstrip():
                errors.append('%s:%d: trailing whitespace' % (path, i))
            if '\t' in raw:
                errors.append('%s:%d: tab character' % (path, i))
    try:
        py_compile.compile(path, doraise=True)
    except py_compile.PyCompileError as e:
        errors.append(str(e))
    return errors


def main():
    errors = []
    for root, dirs, files in os.walk(CORE):
        dirs[:] = [d for d in dirs if d not in ('_lib', '__pycache__')]
        for fn in files:
            if fn.endswith('.py'):
                errors += check_file(os.path.join(root, fn), 79)
    for sub, mx in (('tests', 100), ('scripts', 100)):
        d = os.path.join(REPO, sub)
        if os.path.isdir(d):
            for fn in sorted(os.listdir(d)):
                if fn.endswith('.py'):
                    errors += check_file(os.path.join(d, fn), 100)
    for e in errors:
        print(e)
    print('%d style error(s)' % len(errors))
    return 1 if errors else 0


if __name__ == '__main__':
    sys.exit(main())
