/* Built-in library self-test (replaces reference src/bifrost/testsuite.h:6).
 * Returns the number of failed checks; 0 means the library is healthy. */
#ifndef BF_TESTSUITE_H_INCLUDE_GUARD_
#define BF_TESTSUITE_H_INCLUDE_GUARD_

#ifdef __cplusplus
extern "C" {
#endif

int bfTestSuite(void);

#ifdef __cplusplus
}
#endif

#endif /* BF_TESTSUITE_H_INCLUDE_GUARD_ */
