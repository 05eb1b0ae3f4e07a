/* mname shim: just the Protocol constant tables the reference's dns
 * tests use to build parsed-packet objects by hand. */
'use strict';

module.exports = {
	Protocol: {
		opCodes: { QUERY: 0, IQUERY: 1, STATUS: 2 },
		rCodes: { NOERROR: 0, FORMERR: 1, SERVFAIL: 2,
		    NXDOMAIN: 3, NOTIMP: 4, REFUSED: 5 },
		queryTypes: { A: 1, NS: 2, CNAME: 5, SOA: 6, PTR: 12,
		    MX: 15, TXT: 16, AAAA: 28, SRV: 33, OPT: 41 },
		qClasses: { IN: 1, CS: 2, CH: 3, HS: 4, ANY: 255 }
	}
};
