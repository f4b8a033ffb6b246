import importlib.util


def package_available(name: str) -> bool:
    try:
        return importlib.util.find_spec(name) is not None
    except (ImportError, ValueError, ModuleNotFoundError):
        return False


class RequirementCache:
    def __init__(self, requirement: str = "", module: str = None):
        self.requirement = requirement
        self.module = module or requirement.split(">")[0].split("=")[0].split("<")[0].strip()

    def __bool__(self):
        return package_available(self.module)

    def __str__(self):
        return f"RequirementCache({self.requirement})"

    __repr__ = __str__
