/* shim: this file shadows the kernel-uapi <linux/errno.h> that glibc's
 * <errno.h> includes via <bits/errno.h>, so it must define the errno
 * constants itself (values = asm-generic/errno-base.h ABI). */
#ifndef ROCNR_SHIM_LINUX_ERRNO_H_
#define ROCNR_SHIM_LINUX_ERRNO_H_

#ifndef EPERM
#define EPERM 1
#endif
#ifndef ENOENT
#define ENOENT 2
#endif
#ifndef EIO
#define EIO 5
#endif
#ifndef E2BIG
#define E2BIG 7
#endif
#ifndef EAGAIN
#define EAGAIN 11
#endif
#ifndef ENOMEM
#define ENOMEM 12
#endif
#ifndef EACCES
#define EACCES 13
#endif
#ifndef EFAULT
#define EFAULT 14
#endif
#ifndef EBUSY
#define EBUSY 16
#endif
#ifndef ENODEV
#define ENODEV 19
#endif
#ifndef EINVAL
#define EINVAL 22
#endif
#ifndef ENOTTY
#define ENOTTY 25
#endif
#ifndef ERANGE
#define ERANGE 34
#endif
#ifndef ENOSYS
#define ENOSYS 38
#endif

#include "../rocnr_shim_all.h"

#endif /* ROCNR_SHIM_LINUX_ERRNO_H_ */
