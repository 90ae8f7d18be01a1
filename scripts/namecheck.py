"""Crude undefined-name detector (pyflakes stand-in; none is installed in
this offline image): flags Name loads not bound in any enclosing scope, not
imported and not builtins.  Exists because GPU-only code paths cannot be
executed in CPU CI — a typo there would only surface on the GPU box.
Run: python scripts/namecheck.py <dir>"""
import ast
import builtins
import pathlib
import sys

BUILTINS = set(dir(builtins)) | {"__file__", "__name__", "__doc__",
                                 "__package__", "__spec__", "__loader__",
                                 "__builtins__"}

class Scope:
    def __init__(self, parent=None):
        self.parent = parent
        self.bound = set()
    def bind(self, name): self.bound.add(name)
    def has(self, name):
        s = self
        while s:
            if name in s.bound: return True
            s = s.parent
        return False

def collect_bindings(node, scope):
    # pre-pass: all assignments/defs in this scope body (python hoists names)
    for child in ast.walk(node):
        if isinstance(child, (ast.FunctionDef, ast.AsyncFunctionDef, ast.ClassDef)):
            if child is not node:
                continue
    return scope

class Checker(ast.NodeVisitor):
    def __init__(self, path):
        self.path = path
        self.problems = []
        self.scopes = [Scope()]
    def scope(self): return self.scopes[-1]
    def bind_targets(self, t):
        if isinstance(t, ast.Name): self.scope().bind(t.id)
        elif isinstance(t, (ast.Tuple, ast.List)):
            for e in t.elts: self.bind_targets(e)
        elif isinstance(t, ast.Starred): self.bind_targets(t.value)
    def prebind(self, body):
        for st in body:
            for n in ast.walk(st):
                if isinstance(n, (ast.FunctionDef, ast.AsyncFunctionDef, ast.ClassDef)):
                    self.scope().bind(n.name)
                elif isinstance(n, ast.Assign):
                    for t in n.targets: self.bind_targets(t)
                elif isinstance(n, (ast.AugAssign, ast.AnnAssign)):
                    self.bind_targets(n.target)
                elif isinstance(n, (ast.For, ast.AsyncFor)):
                    self.bind_targets(n.target)
                elif isinstance(n, (ast.Import, ast.ImportFrom)):
                    for a in n.names:
                        self.scope().bind((a.asname or a.name).split('.')[0])
                elif isinstance(n, ast.With):
                    for item in n.items:
                        if item.optional_vars: self.bind_targets(item.optional_vars)
                elif isinstance(n, ast.ExceptHandler):
                    if n.name: self.scope().bind(n.name)
                elif isinstance(n, (ast.comprehension,)):
                    self.bind_targets(n.target)
                elif isinstance(n, ast.NamedExpr):
                    self.bind_targets(n.target)
                elif isinstance(n, (ast.Global, ast.Nonlocal)):
                    for nm in n.names: self.scope().bind(nm)
    def visit_Module(self, node):
        self.prebind(node.body)
        self.generic_visit(node)
    def _visit_func(self, node):
        self.scope().bind(node.name)
        for d in node.decorator_list: self.visit(d)
        sc = Scope(self.scope())
        self.scopes.append(sc)
        a = node.args
        for arg in a.posonlyargs + a.args + a.kwonlyargs:
            sc.bind(arg.arg)
        if a.vararg: sc.bind(a.vararg.arg)
        if a.kwarg: sc.bind(a.kwarg.arg)
        self.prebind(node.body)
        for st in node.body: self.visit(st)
        self.scopes.pop()
    visit_FunctionDef = _visit_func
    visit_AsyncFunctionDef = _visit_func
    def visit_ClassDef(self, node):
        self.scope().bind(node.name)
        for d in node.decorator_list: self.visit(d)
        for b in node.bases: self.visit(b)
        sc = Scope(self.scopes[0])  # class body sees module scope (approx)
        # also allow names from enclosing function scopes (approximation: all)
        sc.parent = self.scope()
        self.scopes.append(sc)
        self.prebind(node.body)
        for st in node.body: self.visit(st)
        self.scopes.pop()
    def visit_Lambda(self, node):
        sc = Scope(self.scope())
        self.scopes.append(sc)
        a = node.args
        for arg in a.posonlyargs + a.args + a.kwonlyargs: sc.bind(arg.arg)
        if a.vararg: sc.bind(a.vararg.arg)
        if a.kwarg: sc.bind(a.kwarg.arg)
        self.visit(node.body)
        self.scopes.pop()
    def _comp(self, node):
        sc = Scope(self.scope())
        self.scopes.append(sc)
        for gen in node.generators:
            self.bind_targets(gen.target)
        self.generic_visit(node)
        self.scopes.pop()
    visit_ListComp = _comp
    visit_SetComp = _comp
    visit_DictComp = _comp
    visit_GeneratorExp = _comp
    def visit_Name(self, node):
        if isinstance(node.ctx, ast.Load):
            if not self.scope().has(node.id) and node.id not in BUILTINS:
                self.problems.append((self.path, node.lineno, node.id))
        else:
            self.scope().bind(node.id)

def check_tree(root):
    problems = []
    for p in sorted(pathlib.Path(root).rglob('*.py')):
        if '__pycache__' in str(p):
            continue
        c = Checker(str(p))
        c.visit(ast.parse(p.read_text()))
        problems.extend(c.problems)
    return problems


if __name__ == '__main__':
    found = check_tree(sys.argv[1])
    for path, line, name in found:
        print(f'{path}:{line}: undefined name {name!r}')
    sys.exit(1 if found else 0)
