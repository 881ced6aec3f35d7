/*
 * Minimal static `rm` for the distroless CC-manager image.
 *
 * The runtime image has no shell and no coreutils, but the image build
 * needs a cleanup step and the DaemonSet's preStop hook must delete the
 * readiness file (same role as the reference's rmsrc/rm.c; this is an
 * independent implementation: recursive descent over openat/fdopendir
 * rather than nftw, so it is immune to path-length limits and does not
 * follow symlinks).
 *
 * Usage: rm [-r] [-f] PATH...
 *   -r  remove directories recursively
 *   -f  ignore missing files and errors
 */

#define _GNU_SOURCE
#include <dirent.h>
#include <errno.h>
#include <fcntl.h>
#include <stdio.h>
#include <string.h>
#include <sys/stat.h>
#include <unistd.h>

static int opt_recursive = 0;
static int opt_force = 0;
static int status = 0;

static void complain(const char *path, const char *what) {
  if (!opt_force) {
    fprintf(stderr, "rm: %s '%s': %s\n", what, path, strerror(errno));
    status = 1;
  }
}

/* Remove everything inside the directory open at dfd, then nothing else
 * (the caller removes the directory entry itself). */
static int empty_dir(int dfd, const char *path) {
  DIR *d = fdopendir(dfd);
  if (!d) {
    close(dfd);
    return -1;
  }
  struct dirent *e;
  int rc = 0;
  while ((e = readdir(d)) != NULL) {
    if (strcmp(e->d_name, ".") == 0 || strcmp(e->d_name, "..") == 0)
      continue;
    if (unlinkat(dirfd(d), e->d_name, 0) == 0)
      continue;
    if (errno != EISDIR && errno != EPERM) {
      complain(e->d_name, "cannot remove");
      rc = -1;
      continue;
    }
    /* descend */
    int sub = openat(dirfd(d), e->d_name, O_RDONLY | O_DIRECTORY | O_NOFOLLOW);
    if (sub < 0) {
      complain(e->d_name, "cannot open");
      rc = -1;
      continue;
    }
    if (empty_dir(sub, e->d_name) != 0)
      rc = -1;
    if (unlinkat(dirfd(d), e->d_name, AT_REMOVEDIR) != 0) {
      complain(e->d_name, "cannot rmdir");
      rc = -1;
    }
  }
  closedir(d); /* also closes dfd */
  (void)path;
  return rc;
}

static void remove_path(const char *path) {
  if (unlink(path) == 0)
    return;
  if (errno == ENOENT) {
    if (!opt_force) {
      fprintf(stderr, "rm: cannot remove '%s': %s\n", path, strerror(errno));
      status = 1;
    }
    return;
  }
  if (errno != EISDIR && errno != EPERM) {
    complain(path, "cannot remove");
    return;
  }
  if (!opt_recursive) {
    errno = EISDIR;
    complain(path, "is a directory");
    return;
  }
  int dfd = open(path, O_RDONLY | O_DIRECTORY | O_NOFOLLOW);
  if (dfd < 0) {
    complain(path, "cannot open");
    return;
  }
  if (empty_dir(dfd, path) != 0 && !opt_force)
    status = 1;
  if (rmdir(path) != 0)
    complain(path, "cannot rmdir");
}

int main(int argc, char **argv) {
  int i = 1;
  for (; i < argc && argv[i][0] == '-' && argv[i][1]; i++) {
    if (strcmp(argv[i], "--") == 0) {
      i++;
      break;
    }
    for (const char *f = argv[i] + 1; *f; f++) {
      switch (*f) {
      case 'r':
      case 'R':
        opt_recursive = 1;
        break;
      case 'f':
        opt_force = 1;
        break;
      default:
        fprintf(stderr, "rm: unknown option -%c\n", *f);
        return 2;
      }
    }
  }
  if (i >= argc) {
    if (opt_force)
      return 0;
    fprintf(stderr, "usage: rm [-rf] PATH...\n");
    return 2;
  }
  for (; i < argc; i++)
    remove_path(argv[i]);
  return status;
}
