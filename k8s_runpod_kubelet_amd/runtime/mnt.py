"""mount(2) flag constants (no python binding exposes them) + helpers for
assembling the child-side mount plan consumed by the native launcher."""

MS_RDONLY = 0x1
MS_NOSUID = 0x2
MS_NODEV = 0x4
MS_NOEXEC = 0x8
MS_BIND = 0x1000
MS_REC = 0x4000
MS_PRIVATE = 0x40000

# (src, dst, fstype, data, flags, readonly) — the native tuple shape
Mount = tuple


def bind(src: str, dst: str, ro: bool = False, rec: bool = True) -> Mount:
    flags = MS_BIND | (MS_REC if rec else 0)
    return (src, dst, "", "", flags, ro)


def proc(dst: str) -> Mount:
    return ("proc", dst, "proc", "", MS_NOSUID | MS_NODEV | MS_NOEXEC, False)


def tmpfs(dst: str, opts: str = "mode=1777") -> Mount:
    return ("tmpfs", dst, "tmpfs", opts, MS_NOSUID | MS_NODEV, False)


def overlay(dst: str, lower: str, upper: str, work: str,
            ro: bool = False) -> Mount:
    data = f"lowerdir={lower},upperdir={upper},workdir={work}"
    return ("overlay", dst, "overlay", data, 0, ro)
